"""Kernel-parameter sweep: run the ablation under each prebuilt variant
(.so dirs under ops/_build_*) in separate processes."""
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.join(HERE, "..")
OPS = os.path.join(ROOT, "dampr_amd", "ops")

variants = [("base", "_build")] + [
    (d[len("_build_"):], d) for d in sorted(os.listdir(OPS))
    if d.startswith("_build_")]

for name, d in variants:
    env = dict(os.environ)
    env["DAMPR_HIP_BUILD_DIR"] = os.path.join(OPS, d)
    print("=== variant", name, flush=True)
    subprocess.run([sys.executable,
                    os.path.join(ROOT, "benchmarks", "ablate_tfidf.py")],
                   env=env, timeout=420)
