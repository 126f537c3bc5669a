"""CPU checks of the C-ABI library: it loads and exports every symbol
include/bydb_gpu.h declares (no compute calls without a GPU)."""
import ctypes
import os
import re

import pytest

import banyandb_amd

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "bydb_gpu.h")
SO = os.path.join(REPO, "banyandb_amd", "libbydb_gpu.so")


def declared_symbols():
    syms = []
    text = open(HEADER).read()
    # function declarations: return type then bydb_... (
    for m in re.finditer(r"\b(bydb_[a-z0-9_]+)\s*\(", text):
        name = m.group(1)
        if name not in syms:
            syms.append(name)
    return syms


def test_so_exports_all_header_symbols():
    lib = ctypes.CDLL(SO)
    missing = []
    for sym in declared_symbols():
        try:
            getattr(lib, sym)
        except AttributeError:
            missing.append(sym)
    assert not missing, f"missing exports: {missing}"


def test_struct_layouts_match_header():
    # ctypes mirrors must track the header structs
    assert ctypes.sizeof(banyandb_amd.BlockDesc) == 144
    assert ctypes.sizeof(banyandb_amd.Partial) == 48
    assert ctypes.sizeof(banyandb_amd.Result) == 72


def test_part_builder_runs_on_cpu():
    b = banyandb_amd.PartBuilder()
    b.gen_series_i64(0, 100, 10 ** 18, 10 ** 6, 0, 1, 7)
    assert b.n_blocks == 1
    assert b.payload_len > 0


def test_session_fails_loudly_without_gpu():
    import torch
    if torch.cuda.is_available():
        return  # covered by GPU tests
    try:
        banyandb_amd.Session(0)
        raise AssertionError("Session() must fail without a GPU")
    except RuntimeError as e:
        assert "no CPU fallback" in str(e) or "failed" in str(e)


def test_header_compiles_as_c99():
    """The drop-in header must stay consumable by cgo: compile and link
    the pure-C consumer (tools/cabi_check.c) with gcc -std=c99."""
    import subprocess, os, tempfile
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    with tempfile.TemporaryDirectory() as td:
        out = os.path.join(td, "cabi_check")
        subprocess.run(
            ["gcc", "-std=c99", "-Wall", "-Werror", "-I",
             os.path.join(repo, "include"),
             os.path.join(repo, "tools", "cabi_check.c"),
             "-L", os.path.join(repo, "banyandb_amd"), "-lbydb_gpu",
             "-Wl,-rpath," + os.path.join(repo, "banyandb_amd"),
             "-o", out],
            check=True)


@pytest.fixture(scope="module")
def cabi_binary(tmp_path_factory):
    import subprocess
    td = tmp_path_factory.mktemp("cabi")
    out = os.path.join(str(td), "cabi_check")
    subprocess.run(
        ["gcc", "-std=c99", "-Wall", "-Werror", "-I",
         os.path.join(REPO, "include"),
         os.path.join(REPO, "tools", "cabi_check.c"),
         "-L", os.path.join(REPO, "banyandb_amd"), "-lbydb_gpu",
         "-Wl,-rpath," + os.path.join(REPO, "banyandb_amd"),
         "-o", out], check=True)
    return out


@pytest.mark.gpu
def test_cabi_consumer_runs_on_gpu(cabi_binary):
    """The pure-C consumer exercises the WHOLE documented surface on a
    real GPU: scalar fold, predicates, per-row group-by, Map partials +
    Combine, frame egress + bydb_reduce_frames dedup, BatchTop, on-disk
    part round trip (the cgo-fidelity check)."""
    import subprocess
    p = subprocess.run([cabi_binary], capture_output=True, text=True,
                       timeout=300)
    print(p.stdout, p.stderr)
    assert p.returncode == 0, p.stderr + p.stdout
    assert "all 7 sections OK" in p.stdout
