#!/bin/bash
# ref_extract.sh — extract the reference's own hot-path scanner code,
# verbatim, from where it lies under /root/reference into
# oracle/_ref/gen/ (gitignored: reference SOURCES never enter this
# repo's history; the committed artifact is this recipe).
#
# Extracted regions (validated by signature greps below):
#   gamma_index_ivfpq.h:76-389   QueryTables (table builders for
#                                use_precomputed_table 0/1/2, L2 + IP)
#                                + WrappedSearchResult
#   gamma_index_ivfpq.h:409-421  IVFPQScannerT::dis0 + init_list
#   gamma_index_ivfpq.h:923-953  the Gamma ADC scan loop
#                                (scan_list_with_table: delete mask,
#                                IsValid, m-ordered table adds)
#   gamma_index_flat.cc:48-130   FlatScanCtx + ComputeScoreBatch +
#                                FlatScanRange (FLAT scoring + heap)
#   gamma_index_ivfflat.h:36-92  GammaIVFFlatScanner (IVFFLAT scan)
set -euo pipefail

REF=${REF:-/root/reference}
OUT=$(dirname "$0")/_ref/gen
IMPL=$REF/internal/engine/index/impl

if [ ! -d "$IMPL" ]; then
  echo "reference not present at $REF — skipping extraction" >&2
  exit 2
fi

mkdir -p "$OUT"

need() { # need <file> <pattern> — refuse to extract from a drifted ref
  grep -q "$2" "$1" || { echo "MISMATCH: $2 not in $1" >&2; exit 1; }
}

H=$IMPL/gamma_index_ivfpq.h
need "$H" "^struct QueryTables {"
need "$H" "void scan_list_with_table(size_t ncode, const uint8_t \*codes,"
sed -n '76,389p'  "$H" > "$OUT/qtables.inc"
sed -n '409,421p' "$H" > "$OUT/ivfpq_init_list.inc"
sed -n '923,953p' "$H" > "$OUT/gamma_scan.inc"
grep -q "struct QueryTables"          "$OUT/qtables.inc"
grep -q "struct WrappedSearchResult"  "$OUT/qtables.inc"
grep -q "precompute_list_tables_L2"   "$OUT/qtables.inc"
grep -q "void init_list"              "$OUT/ivfpq_init_list.inc"
grep -q "scan_list_with_table"        "$OUT/gamma_scan.inc"
grep -q "kDelIdxMask"                 "$OUT/gamma_scan.inc"
# the extracted regions must be brace-balanced translation units
for f in qtables.inc gamma_scan.inc; do
  python3 - "$OUT/$f" <<'EOF'
import sys
s = open(sys.argv[1]).read()
assert s.count('{') == s.count('}'), f"unbalanced braces in {sys.argv[1]}"
EOF
done

F=$IMPL/gamma_index_flat.cc
need "$F" "^struct FlatScanCtx {"
need "$F" "inline void ComputeScoreBatch"
sed -n '48,130p' "$F" > "$OUT/flat_scan.inc"
grep -q "struct FlatScanCtx"   "$OUT/flat_scan.inc"
grep -q "ComputeScoreBatch"    "$OUT/flat_scan.inc"
grep -q "FlatScanRange"        "$OUT/flat_scan.inc"

V=$IMPL/gamma_index_ivfflat.h
need "$V" "struct GammaIVFFlatScanner : faiss::InvertedListScanner {"
sed -n '35,92p' "$V" > "$OUT/ivfflat_scanner.inc"
grep -q "^template" "$OUT/ivfflat_scanner.inc"
grep -q "GammaIVFFlatScanner"  "$OUT/ivfflat_scanner.inc"
grep -q "scan_codes"           "$OUT/ivfflat_scanner.inc"

echo "extracted $(wc -l "$OUT"/*.inc | tail -1 | awk '{print $1}') reference lines into $OUT" >&2
