"""API-compatibility oracle for HorovodRunner.

Asserts the exact contracts the reference test suite locks
(reference tests/horovod/runner_base_test.py:24-59): byte-for-byte
FullArgSpec of __init__/run, keyword-only enforcement, in-process
execution for np=-1, and return-value passthrough.
"""

from inspect import getfullargspec, FullArgSpec
import unittest

from sparkdl import HorovodRunner


class HorovodRunnerBaseTestCase(unittest.TestCase):

    def test_func_signature(self):
        """__init__ and run signatures match the reference oracle."""
        init_spec = getfullargspec(HorovodRunner.__init__)
        self.assertEqual(init_spec, FullArgSpec(
            args=['self'], varargs=None, varkw=None, defaults=None,
            kwonlyargs=['np', 'driver_log_verbosity'],
            kwonlydefaults={'driver_log_verbosity': 'log_callback_only'},
            annotations={}))
        run_spec = getfullargspec(HorovodRunner.run)
        self.assertEqual(run_spec, FullArgSpec(
            args=['self', 'main'], varargs=None, varkw='kwargs',
            defaults=None, kwonlyargs=[], kwonlydefaults=None,
            annotations={}))

    def test_init_keyword_only(self):
        """np must be passed as a keyword argument."""
        with self.assertRaises(TypeError):
            HorovodRunner(2)

    def test_run(self):
        """np=-1 invokes main in the same process (side effect visible)."""
        hr = HorovodRunner(np=-1)
        data = []

        def append(value):
            data.append(value)

        hr.run(append, value=1)
        self.assertEqual(data[0], 1)

    def test_return_value(self):
        """The return value is returned to the user."""
        hr = HorovodRunner(np=-1)
        return_value = hr.run(lambda: 42)
        self.assertEqual(return_value, 42)

    def test_version(self):
        import sparkdl
        self.assertEqual(sparkdl.__version__, '2.2.0-db1')

    def test_bad_verbosity(self):
        with self.assertRaises(ValueError):
            HorovodRunner(np=-1, driver_log_verbosity="everything")


if __name__ == "__main__":
    unittest.main()
