"""Build: `python setup.py build_ext --inplace` compiles the gfx950 HIP
extension in-tree (equivalent to __graft_entry__.build())."""
import os

from setuptools import find_packages, setup


class _BuildExtCommand:
    pass


def _build_inplace():
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from sparkflow_amd.ops import loader

    loader.build(verbose=True)


if __name__ == "__main__":
    import sys

    if "build_ext" in sys.argv:
        _build_inplace()
        sys.argv = [a for a in sys.argv if a not in ("build_ext", "--inplace")]
        if len(sys.argv) == 1:
            sys.exit(0)

    setup(
        name="sparkflow-amd",
        version="0.1.0",
        description=("MI355X-native asynchronous parameter-server training "
                     "framework with SparkFlow's Spark-ML API"),
        packages=find_packages(include=["sparkflow_amd", "sparkflow_amd.*"]),
        python_requires=">=3.8",
        install_requires=["numpy", "dill"],
        extras_require={"spark": ["pyspark>=2.4"]},
    )
