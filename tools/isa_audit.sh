#!/bin/bash
# Static ISA audit of a HIP kernel (no GPU needed): compiles the source
# to gfx950 assembly and histograms instruction classes for a named
# kernel symbol.  Used to diagnose issue-bound kernels (see
# profiles/r01_chunk_kernel_notes.md §Static ISA audit).
#
# usage: tools/isa_audit.sh <file.hip> <symbol-substring>
set -e
SRC=${1:?usage: isa_audit.sh <file.hip> <symbol-substring>}
SYM=${2:?kernel symbol substring}
HIPDIR=$(dirname "$SRC")
OUT=/tmp/isa_audit_$$.s
TORCH_INC=$(python3 -c "import torch.utils.cpp_extension as c; print(' '.join('-I'+p for p in c.include_paths()))")
/opt/rocm/bin/hipcc --offload-arch=gfx950 -O3 -S --offload-device-only \
  "$SRC" -o "$OUT" -I"$HIPDIR" -I/usr/include/python3.10 $TORCH_INC \
  -D__HIP_PLATFORM_AMD__ 2>/dev/null
LABEL=$(grep -oE "^_ZN[0-9a-zA-Z_]*${SYM}[0-9a-zA-Z_]*:" "$OUT" | head -1 | tr -d ':')
if [ -z "$LABEL" ]; then echo "symbol not found: $SYM"; exit 1; fi
START=$(grep -n "^${LABEL}:" "$OUT" | head -1 | cut -d: -f1)
BODY=$(awk -v s="$START" 'NR>=s && /s_endpgm/{exit} NR>=s' "$OUT")
TOTAL=$(echo "$BODY" | grep -cE "^\s+[a-z]")
echo "kernel: $LABEL"
echo "static instructions: $TOTAL"
echo "$BODY" | grep -oE "^\s+[a-z][a-z0-9_]+" | sort | uniq -c | sort -rn | head -25
rm -f "$OUT"
