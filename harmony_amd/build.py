"""Build libhbls.so (in-tree, so the .so travels with the repo snapshot).

  python -m harmony_amd.build

hipcc cross-compiles for gfx950 without a GPU present."""
import os
import subprocess
import sys

DIR = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(DIR, "csrc", "hbls_dev.hip")
OUT = os.path.join(DIR, "libhbls.so")

CMD = [
    "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-w",
    "-shared", "-fPIC", SRC, "-o", OUT,
]
# instrumented twin (fp_mul call counter; loaded explicitly by experiments)
OUT_COUNT = os.path.join(DIR, "libhbls_count.so")
CMD_COUNT = CMD[:-1] + [OUT_COUNT, "-DHBLS_COUNT_MULS"]
# compact-code twin: fp2 ops call the register-ABI CIOS instead of inlining it
OUT_CS = os.path.join(DIR, "libhbls_cs.so")
CMD_CS = CMD[:-1] + [OUT_CS, "-DHBLS_FP2_RS"]


def build_cs(force=False):
    if not force and os.path.exists(OUT_CS) and \
            os.path.getmtime(OUT_CS) > os.path.getmtime(SRC):
        return OUT_CS
    print("+", " ".join(CMD_CS), flush=True)
    subprocess.check_call(CMD_CS)
    return OUT_CS


def build_count(force=False):
    if not force and os.path.exists(OUT_COUNT) and \
            os.path.getmtime(OUT_COUNT) > os.path.getmtime(SRC):
        return OUT_COUNT
    print("+", " ".join(CMD_COUNT), flush=True)
    subprocess.check_call(CMD_COUNT)
    return OUT_COUNT


def build(force=False):
    if not force and os.path.exists(OUT) and \
            os.path.getmtime(OUT) > os.path.getmtime(SRC):
        return OUT
    print("+", " ".join(CMD), flush=True)
    subprocess.check_call(CMD)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print("built", OUT)
