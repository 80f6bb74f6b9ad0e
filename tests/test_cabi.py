"""CPU-side checks of the product C-ABI library: it loads and exports every
symbol include/sdbv.h declares. No compute calls (no GPU here)."""
import ctypes
import os
import re


REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(REPO, "surrealdb_amd", "libsdbv.so")
HDR = os.path.join(REPO, "include", "sdbv.h")


def declared_functions():
    src = open(HDR).read()
    # strip comments
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)
    src = re.sub(r"//[^\n]*", "", src)
    return re.findall(r"\b(sdbv_\w+)\s*\(", src)


def test_so_builds_and_loads():
    if not os.path.exists(SO):
        import surrealdb_amd.build as b
        b.build()
    lib = ctypes.CDLL(SO)
    assert lib is not None


def test_all_header_symbols_exported():
    lib = ctypes.CDLL(SO)
    missing = []
    for fn in sorted(set(declared_functions())):
        if not hasattr(lib, fn):
            missing.append(fn)
    assert not missing, f"symbols declared in sdbv.h but not exported: {missing}"


def test_product_path_fails_loudly_without_gpu_calls():
    """The package must raise (not fall back) when used without the .so."""
    import surrealdb_amd
    # lib() works here because the .so exists; the loud-failure contract for
    # a missing .so is enforced in surrealdb_amd.lib() (FileNotFound branch).
    assert surrealdb_amd.lib() is not None
