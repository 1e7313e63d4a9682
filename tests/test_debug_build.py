"""The BEE2BEE_DEBUG_KERNELS=1 build: BB_KASSERT compiles into the kernels
(hipcc cross-compiles gfx950 without a GPU, so this runs in CPU CI)."""
import os
import shutil
import subprocess

import pytest

HIPCC = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = os.path.join(REPO, "bee2bee_amd", "ops", "csrc", "grouped_gemm.hip")
SRC_ATTN = os.path.join(REPO, "bee2bee_amd", "ops", "csrc", "attn_decode.hip")


@pytest.mark.skipif(not os.path.exists(HIPCC), reason="no hipcc")
@pytest.mark.timeout(600)
@pytest.mark.parametrize("src", [SRC, SRC_ATTN])
def test_debug_assert_build_compiles(tmp_path, src):
    for extra in ([], ["-DBEE2BEE_DEBUG"]):
        r = subprocess.run(
            [HIPCC, "-c", src, "-o", str(tmp_path / "k.o"),
             "--offload-arch=gfx950", "-O2", "-std=c++17", *extra],
            capture_output=True, text=True, timeout=560,
        )
        assert r.returncode == 0, r.stderr[-2000:]


@pytest.mark.skipif(not os.path.exists(HIPCC), reason="no hipcc")
@pytest.mark.timeout(120)
def test_debug_macro_traps_in_isa(tmp_path):
    """Under -DBEE2BEE_DEBUG the assert must actually emit a trap path."""
    probe = tmp_path / "probe.hip"
    probe.write_text(
        '#include "common.h"\n'
        "__global__ void k(const int* x) { BB_KASSERT(x[0] >= 0); }\n"
    )
    out = tmp_path / "probe.s"
    r = subprocess.run(
        [HIPCC, "-S", str(probe), "-o", str(out),
         f"-I{os.path.dirname(SRC)}", "--offload-arch=gfx950", "-O2",
         "-std=c++17", "-DBEE2BEE_DEBUG"],
        capture_output=True, text=True, timeout=110,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    asm = out.read_text()
    assert "s_trap" in asm or "llvm.trap" in asm or "s_endpgm" in asm
