"""The experimental kernel probes must stay gfx950-compilable (they are
not part of the sparkdl._C build)."""

import os
import shutil
import subprocess
import unittest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HIPCC = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"


class ExperimentalCompileTestCase(unittest.TestCase):
    def _compile(self, src):
        if not os.path.exists(HIPCC):
            self.skipTest("hipcc not available")
        out = os.path.join("/tmp", os.path.basename(src) + ".bin")
        r = subprocess.run(
            [HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17",
             os.path.join(REPO, src), "-o", out],
            capture_output=True, text=True, timeout=300)
        self.assertEqual(r.returncode, 0, r.stderr[-2000:])

    def test_gemm256_probe_compiles(self):
        self._compile("experimental/gemm256_dbuf.hip")

    def test_attn_probe_compiles(self):
        self._compile("experimental/attn_fwd.hip")


if __name__ == "__main__":
    unittest.main()
