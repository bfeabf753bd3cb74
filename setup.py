"""Install dampr_amd (API parity with the reference's setup.py; the
reference installs a pure-Python package, this one additionally carries
gfx950 HIP sources).

The HIP extension builds IN-TREE on first use (dampr_amd/ops/native.py)
or explicitly via:

    python -c "import __graft_entry__ as g; g.build()"

so the compiled .so lives under dampr_amd/ops/_build and travels with the
source tree; setup.py deliberately does not relocate it into
site-packages.
"""
from setuptools import find_packages, setup

setup(
    name="dampr_amd",
    version="0.1.0",
    description="MI355X-native out-of-core dataflow engine with the "
                "Dampr API",
    packages=find_packages(include=["dampr_amd", "dampr_amd.*"]),
    package_data={"dampr_amd.ops": ["hip/*.hip", "hip/*.h"]},
    python_requires=">=3.8",
    test_suite="tests",
)
