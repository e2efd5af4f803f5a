#!/usr/bin/env bash
# Cross-check tests/golden/oracle_kats.json against the REAL harmony-one
# libbls (herumi bls + mcl forks, cgo), the library the reference node links
# (reference Makefile:71-73 `make -C bls BLS_SWAP_G=1`, Dockerfile:34-40).
#
# This is the one-command closure of the round-1 parity gap (SURVEY.md §8c,
# DESIGN.md §2): the reference ships no sig/hash KATs, so the Fp2 sqrt root
# choice and the default G2 cofactor-clearing mode of the harmony-one/mcl
# fork are pinned only by restatement.  Run this the moment a network + Go
# toolchain exist; it reports, per KAT category and per cofactor mode,
# whether the committed vectors are bit-exact against the real cgo path.
#
# Requirements (NOT present in the build container — that is the point):
#   go >= 1.16, gcc/g++, make, git, network access (or pre-cloned repos).
#
# Usage:
#   tools/crosscheck_libbls.sh [WORKDIR]
# Environment:
#   BLS_REPO / MCL_REPO   override clone URLs (or point at local clones)
#   BLS_REF               git ref for harmony-one/bls (default v0.0.6, the
#                         go.mod pin — reference go.mod:27)
#   MCL_REF               git ref for harmony-one/mcl (default master: the
#                         reference Dockerfile clones master, unpinned)
set -euo pipefail

HERE="$(cd "$(dirname "$0")" && pwd)"
REPO_ROOT="$(dirname "$HERE")"
KATS="$REPO_ROOT/tests/golden/oracle_kats.json"
GOLDEN_SK_PK="$REPO_ROOT/tests/golden/sk_pk.json"
WORK="${1:-$REPO_ROOT/gpurun_out/crosscheck}"
BLS_REPO="${BLS_REPO:-https://github.com/harmony-one/bls.git}"
MCL_REPO="${MCL_REPO:-https://github.com/harmony-one/mcl.git}"
BLS_REF="${BLS_REF:-v0.0.6}"
MCL_REF="${MCL_REF:-master}"

[ -f "$KATS" ] || { echo "missing $KATS" >&2; exit 1; }
command -v go >/dev/null || { echo "go toolchain required" >&2; exit 1; }

mkdir -p "$WORK"
cd "$WORK"

# 1. clone + pin
[ -d mcl ] || git clone "$MCL_REPO" mcl
[ -d bls ] || git clone "$BLS_REPO" bls
git -C mcl checkout -q "$MCL_REF"
git -C bls checkout -q "$BLS_REF"

# 2. build exactly like the reference (Makefile `libs` target)
make -C mcl -j"$(nproc)"
make -C bls BLS_SWAP_G=1 -j"$(nproc)"

# 3. build + run the Go diff program against the committed KATs
mkdir -p gocheck && cp "$HERE/crosscheck_libbls.go" gocheck/main.go
cd gocheck
[ -f go.mod ] || {
    go mod init crosscheck
    go mod edit -require=github.com/harmony-one/bls@"$BLS_REF"
    go mod edit -replace=github.com/harmony-one/bls="$WORK/bls"
    go mod tidy
}
export CGO_CFLAGS="-I$WORK/bls/include -I$WORK/mcl/include"
export CGO_LDFLAGS="-L$WORK/bls/lib -L$WORK/mcl/lib"
export LD_LIBRARY_PATH="$WORK/bls/lib:$WORK/mcl/lib:${LD_LIBRARY_PATH:-}"
go build -o crosscheck .
./crosscheck -kats "$KATS" ${GOLDEN_SK_PK:+-skpk "$GOLDEN_SK_PK"} | tee "$WORK/crosscheck_report.txt"

# Expected-diff format (what the report prints per category):
#   sk_pk:       OK n=…            every committed sk->pk matches the cgo path
#   hash_to_g2:  OK mode=fast|full  which committed cofactor-mode column the
#                                   real fork's SignHash(sk=1) output equals —
#                                   this single line settles the G2 cofactor
#                                   default AND the Fp2 sqrt convention
#   signatures:  OK n=…            SignHash bit-exact + VerifyHash accepts
#   aggregate:   OK                aggregate sig + masked-key verify agrees
# Any MISMATCH line prints the index, the committed hex and the cgo hex.
# If hash_to_g2 matches the `full_h2` column instead of `fast`, flip the
# default with hbls_set_g2_cofactor_mode(0) / oracle_set_g2_cofactor_mode(0)
# and re-run the repo's parity suite: both modes are implemented and tested.
