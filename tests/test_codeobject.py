"""Static checks on the SHIPPED gfx950 code object (CPU-side, no GPU).

Extracts the gfx950 hsaco from the in-tree extension's fat binary and
disassembles it: the GEMM kernels must actually be MFMA kernels (matrix
instructions present, LDS-staged, barriered) — a guard against any silent
fallback to scalar code, complementing the GPU-side numerics tests.
"""

import os
import re
import struct
import subprocess

import pytest

SO_PATH = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                       "sparse_coding_amd", "ops", "_sae_hip.so")
OBJDUMP = "/opt/rocm/lib/llvm/bin/llvm-objdump"


def _extract_hsaco(tmp_path):
    data = open(SO_PATH, "rb").read()
    magic = b"__CLANG_OFFLOAD_BUNDLE__"
    i = data.find(magic)
    assert i >= 0, "no offload bundle in _sae_hip.so"
    off = i + len(magic)
    (num,) = struct.unpack_from("<Q", data, off)
    off += 8
    for _ in range(num):
        o, sz, idl = struct.unpack_from("<QQQ", data, off)
        off += 24
        ident = data[off:off + idl].decode()
        off += idl
        if "gfx950" in ident and sz > 0:
            p = tmp_path / "k950.hsaco"
            p.write_bytes(data[i + o:i + o + sz])
            return str(p)
    raise AssertionError("no gfx950 code object found in the fat binary")


@pytest.fixture(scope="module")
def disasm(tmp_path_factory):
    if not os.path.exists(SO_PATH):
        pytest.skip("extension not built (run python -m sparse_coding_amd.ops.build)")
    if not os.path.exists(OBJDUMP):
        pytest.skip("llvm-objdump not available")
    hsaco = _extract_hsaco(tmp_path_factory.mktemp("cobj"))
    out = subprocess.run([OBJDUMP, "-d", "--mcpu=gfx950", hsaco],
                         capture_output=True, text=True, check=True).stdout
    # split per kernel symbol
    sections = {}
    cur = None
    for line in out.splitlines():
        m = re.match(r"^[0-9a-f]+ <(.+)>:$", line)
        if m:
            cur = m.group(1)
            sections[cur] = []
        elif cur is not None:
            sections[cur].append(line)
    return {k: "\n".join(v) for k, v in sections.items()}


GEMM_KERNELS = ["k_enc_fwd_t", "k_dec_fwd_t", "k_gc_t", "k_grad_w_t",
                "k_enc_fwd2_t", "k_gc2_t", "k_gc_thresh_t"]


def _find(disasm, stem):
    return [body for name, body in disasm.items() if stem in name]


@pytest.mark.parametrize("stem", GEMM_KERNELS)
def test_gemm_kernels_are_mfma(disasm, stem):
    bodies = _find(disasm, stem)
    assert len(bodies) >= 2, f"{stem}: both TBK variants must be compiled"
    for body in bodies:
        assert "v_mfma_f32_32x32x2" in body, f"{stem}: no f32 MFMA instructions"
        assert "ds_write" in body and "ds_read" in body, f"{stem}: no LDS staging"
        assert "s_barrier" in body, f"{stem}: no barrier (pipelined staging gone?)"


def test_mfma_count_matches_tiling(disasm):
    """Each K-step of the 128x128 tile issues TBK/2 * 2 MFMA per wave; the
    unrolled loop body must contain at least TBK MFMA instructions."""
    for name, body in disasm.items():
        if "k_enc_fwd_tILi16" in name:
            assert body.count("v_mfma_f32_32x32x2") >= 16
        if "k_enc_fwd_tILi32" in name:
            assert body.count("v_mfma_f32_32x32x2") >= 32


def test_small_kernels_present(disasm):
    for stem in ("k_row_norms", "k_project_adam", "k_bias_adam",
                 "k_transpose_scale", "k_lista_bwd_elem", "k_resample",
                 "k_topk_select"):
        assert _find(disasm, stem), f"{stem} missing from code object"
