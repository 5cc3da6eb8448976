"""Smoke tests: the CPU-runnable examples execute end-to-end."""

import os
import subprocess
import sys
import unittest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class ExamplesTestCase(unittest.TestCase):
    def _run(self, *argv):
        return subprocess.run(
            [sys.executable] + list(argv), cwd=REPO, timeout=600,
            capture_output=True, text=True)

    def test_mnist_mlp_example(self):
        r = self._run("examples/mnist_mlp.py", "--np", "-2",
                      "--epochs", "1")
        self.assertEqual(r.returncode, 0, r.stdout + r.stderr)
        self.assertIn("rank-0 returned final loss", r.stdout)
        self.assertIn("Epoch 0", r.stdout)  # LogCallback via log_to_driver

    def test_xgboost_example(self):
        r = self._run("examples/xgboost_tabular.py")
        self.assertEqual(r.returncode, 0, r.stdout + r.stderr)
        self.assertIn("save/load roundtrip ok", r.stdout)


if __name__ == "__main__":
    unittest.main()
