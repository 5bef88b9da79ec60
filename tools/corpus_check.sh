#!/bin/bash
# corpus_check.sh — one-command external parity pin (INTEGRATION.md §
# "Verifying against real Ceph").
#
# Usage:
#   tools/corpus_check.sh <corpus-dir> [plugin]
#
# <corpus-dir> holds directories in ceph-erasure-code-corpus layout
# (`plugin=X stripe-width=W k=K m=M ...` with chunk files 0..n-1 and
# `content`), e.g. a checkout of ceph/ceph-erasure-code-corpus or a
# directory produced on a real Ceph install with
# ceph_erasure_code_non_regression --create. Every directory whose
# parameters this backend supports is replayed against [plugin]
# (default mi355x — needs a GPU; use `oracle` for the CPU restatement).
#
# Exit 0 = every supported directory re-encodes and re-decodes
# byte-identically; any mismatch prints the failing directory and chunk.
set -u
ROOT="$(cd "$(dirname "$0")/.." && pwd)"
TOOL="$ROOT/ceph_amd/harness/ec_non_regression"
PLUGDIR="$ROOT/ceph_amd/harness"
BASE="${1:?usage: corpus_check.sh <corpus-dir> [plugin]}"
PLUGIN="${2:-mi355x}"

[ -x "$TOOL" ] || { echo "build first: make harness"; exit 2; }

# plugin-name mapping: upstream corpora are written by jerasure/isa/lrc/
# shec/clay plugins; this backend serves jerasure+isa techniques through
# the single 'mi355x' plugin (oracle for CPU checks), and lrc/shec/clay
# through its composite plugins.
map_plugin() {
  case "$1" in
    jerasure|isa) echo "$PLUGIN" ;;
    lrc|shec|clay) echo "$1" ;;
    mi355x|oracle) echo "$PLUGIN" ;;
    *) echo "" ;;
  esac
}

pass=0; fail=0; skip=0
shopt -s nullglob
for dir in "$BASE"/plugin=*; do
  name="$(basename "$dir")"
  src_plugin="${name#plugin=}"; src_plugin="${src_plugin%% *}"
  dst_plugin="$(map_plugin "$src_plugin")"
  if [ -z "$dst_plugin" ]; then
    echo "SKIP $name (no equivalent plugin)"; skip=$((skip+1)); continue
  fi
  args=()
  for tok in $name; do
    case "$tok" in
      plugin=*) ;;
      stripe-width=*) args+=(-s "${tok#stripe-width=}") ;;
      technique=reed_sol_van)
        # name collision: plugin=jerasure's reed_sol_van is the jerasure
        # Vandermonde matrix, served here as jerasure_reed_sol_van; the
        # isa plugin's reed_sol_van keeps its name
        if [ "$src_plugin" = jerasure ]; then
          args+=(-P technique=jerasure_reed_sol_van)
        else
          args+=(-P "$tok")
        fi ;;
      *) args+=(-P "$tok") ;;
    esac
  done
  # the tool keys the directory name on the writing plugin: stage a
  # renamed copy so mi355x reads corpora written by jerasure/isa
  workdir="$BASE"
  if [ "$src_plugin" != "$dst_plugin" ]; then
    workdir="$(mktemp -d)"
    cp -r "$dir" "$workdir/$(echo "$name" | sed "s/^plugin=$src_plugin/plugin=$dst_plugin/")"
  fi
  if "$TOOL" -d "$PLUGDIR" --base "$workdir" -p "$dst_plugin" "${args[@]}" --check; then
    echo "OK   $name"
    pass=$((pass+1))
  else
    echo "FAIL $name"
    fail=$((fail+1))
  fi
  [ "$workdir" != "$BASE" ] && rm -rf "$workdir"
done
echo "corpus check: $pass ok, $fail failed, $skip skipped"
[ "$fail" -eq 0 ]
