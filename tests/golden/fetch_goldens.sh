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
chmod -R u+w "$DST"/legacy_oa_*
