#!/usr/bin/env bash
# Stage mlsl_amd into an install prefix (reference scripts/install.sh
# analog): lib/, include/mlsl/, python package, samples and env script.
set -euo pipefail
PREFIX="${1:-/opt/mlsl_amd}"
ROOT="$(cd "$(dirname "$0")/.." && pwd)"

make -C "$ROOT" -j lib samples apitest

mkdir -p "$PREFIX"/{lib,include/mlsl,python/mlsl_amd,bin,doc}
cp "$ROOT"/mlsl_amd/libmlsl_amd.so            "$PREFIX"/lib/
cp "$ROOT"/mlsl_amd/csrc/include/mlsl/*.h*    "$PREFIX"/include/mlsl/
cp -r "$ROOT"/mlsl_amd/*.py "$ROOT"/mlsl_amd/ops "$ROOT"/mlsl_amd/parallel \
      "$ROOT"/mlsl_amd/models "$ROOT"/mlsl_amd/utils "$PREFIX"/python/mlsl_amd/
cp "$PREFIX"/lib/libmlsl_amd.so               "$PREFIX"/python/mlsl_amd/
cp "$ROOT"/build/mlsl_sample "$ROOT"/build/cmlsl_sample "$ROOT"/build/api_selftest \
      "$PREFIX"/bin/ 2>/dev/null || true
cp "$ROOT"/docs/*.md "$PREFIX"/doc/
sed "s|@PREFIX@|$PREFIX|g" "$ROOT"/scripts/mlslvars.sh > "$PREFIX"/mlslvars.sh

echo "installed to $PREFIX — source $PREFIX/mlslvars.sh to use"
