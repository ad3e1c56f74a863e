"""Invoke-style task entry points (parity with the reference's tasks.py). Usable
without the `invoke` package: `python tasks.py <test|gpu-test|build|bench|check>`."""
import subprocess
import sys


def test():
    return subprocess.call([sys.executable, "-m", "pytest", "tests/", "-x", "-q", "-m", "not gpu"])


def gpu_test():
    return subprocess.call([sys.executable, "-m", "pytest", "tests/", "-x", "-q", "-m", "gpu"])


def build():
    return subprocess.call([sys.executable, "-m", "perceiver_amd.ops.build"])


def bench():
    return subprocess.call([sys.executable, "bench.py", "--steps", "10", "--warmup", "3"])


def check():
    return subprocess.call([sys.executable, "-m", "compileall", "-q", "perceiver_amd", "tests"])


TASKS = {"test": test, "gpu-test": gpu_test, "build": build, "bench": bench, "check": check}

if __name__ == "__main__":
    name = sys.argv[1] if len(sys.argv) > 1 else "test"
    sys.exit(TASKS[name]())
