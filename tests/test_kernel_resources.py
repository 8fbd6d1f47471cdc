"""Static resource audit of the built HIP extension: no kernel may spill to
scratch.

Register spills cost this project twice before they were caught: the
"register-cached" scan kernels spilled at behaviour-batch row counts
(94-148 us on [16384, 1024] LayerNorm backward), and the vectorized and
channels-last kernels spilled 68-342 scratch ops because without
``__launch_bounds__`` the compiler budgets VGPRs for 1024-thread blocks
(128/wave).  llvm-objdump on the gfx950 code object catches both classes
without a GPU, so this runs in the CPU suite.
"""

import re
import shutil
import subprocess
from pathlib import Path

import pytest

LLVM = "/opt/rocm/lib/llvm/bin"
TARGET = "hipv4-amdgcn-amd-amdhsa--gfx950"


def _find_ext():
    ops_dir = Path(__file__).resolve().parent.parent / "sheeprl_amd" / "ops"
    sos = list(ops_dir.glob("_sheep_hip*.so"))
    return sos[0] if sos else None


def test_no_kernel_spills_to_scratch(tmp_path):
    so = _find_ext()
    if so is None:
        pytest.skip("extension not built in-tree")
    if not (Path(LLVM) / "clang-offload-bundler").exists():
        pytest.skip("ROCm LLVM tools unavailable")
    data = so.read_bytes()
    idx = data.find(b"__CLANG_OFFLOAD_BUNDLE__")
    assert idx >= 0, "no offload bundle in the extension"
    bundle = tmp_path / "bundle.bin"
    bundle.write_bytes(data[idx:])
    hsaco = tmp_path / "gfx950.hsaco"
    subprocess.run(
        [f"{LLVM}/clang-offload-bundler", "--unbundle", "--type=o",
         f"--input={bundle}", f"--targets={TARGET}", f"--output={hsaco}"],
        check=True,
    )
    dis = subprocess.run(
        [f"{LLVM}/llvm-objdump", "-d", "--mcpu=gfx950", str(hsaco)],
        check=True, capture_output=True, text=True,
    ).stdout
    cur, count, spilling = None, 0, {}
    for ln in dis.splitlines():
        m = re.match(r"^[0-9a-f]+ <(.+)>:", ln)
        if m:
            if cur and count:
                spilling[cur] = count
            cur, count = m.group(1), 0
        elif "scratch_" in ln:
            count += 1
    if cur and count:
        spilling[cur] = count
    assert not spilling, (
        f"{len(spilling)} kernels use scratch (register spill): "
        + "; ".join(f"{n[:80]}={c}" for n, c in list(spilling.items())[:5])
    )
