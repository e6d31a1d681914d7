#!/bin/bash
# Copies the reference's committed oa-format sstable fixtures into tests/golden/
# so parity tests can run where /root/reference is absent (the GPU box).
# Provenance: apache/cassandra test/data/legacy-sstables/oa/legacy_tables/*,
# committed binary test DATA (not source code), read-verified by the reference's
# own io/sstable/LegacySSTableTest.java.
set -e
SRC=/root/reference/test/data/legacy-sstables/oa/legacy_tables
DST=$(dirname "$0")
for t in legacy_oa_simple legacy_oa_clust; do
  mkdir -p "$DST/$t"
  cp "$SRC/$t"/oa-1-big-* "$DST/$t/"
done
# BTI (version da) fixtures, same provenance (read-verified by the
# reference's LegacySSTableTest): anchors for the round-2 BTI reader/writer
SRCD=/root/reference/test/data/legacy-sstables/da/legacy_tables
for t in legacy_da_simple legacy_da_clust; do
  mkdir -p "$DST/$t"
  cp "$SRCD/$t"/da-1-bti-* "$DST/$t/"
done
chmod -R u+w "$DST"/legacy_oa_* "$DST"/legacy_da_*
# counter-table fixtures (same provenance): anchors for round-2 counter support
for t in legacy_oa_simple_counter legacy_oa_clust_counter; do
  mkdir -p "$DST/$t"
  cp "$SRC/$t"/oa-1-big-* "$DST/$t/"
done
