"""Manual GPU check (round-2 queue): GPU snappy generate parity vs the
oracle generator, then compact+verify of the result. Not yet in the pytest
suite: the last GPU minutes of round 1 ran out before it could be validated
(the writer path it exercises is the same snappy_out writer the green
test_snappy_pipeline already covers)."""
import os, subprocess, sys, tempfile
sys.path.insert(0, os.getcwd())
import cassandra_amd as ca
O = os.path.join(os.getcwd(), "oracle", "bin", "oracle_tool")
CS = ["Data.db","Index.db","CompressionInfo.db","Filter.db","Digest.crc32","Statistics.db","Summary.db","TOC.txt"]
d = tempfile.mkdtemp()
os.makedirs(d+"/o"); os.makedirs(d+"/g")
subprocess.run([O,"gen",d+"/o","seed=11","n=2","rows=2000","vlen=300","overlap=15","tomb=10","snappy=1"],check=True,capture_output=True)
ca.generate(d+"/g", seed=11, n_sstables=2, rows_per_sstable=2000, value_len=300, overlap_pct=15, tombstone_pct=10, snappy=True)
for g in (1,2):
    for c in CS:
        a=open(f"{d}/o/oa-{g}-big-{c}","rb").read(); b=open(f"{d}/g/oa-{g}-big-{c}","rb").read()
        assert a==b, f"gen mismatch oa-{g} {c}"
out=f"{d}/g/oa-90-big"
ca.compact([f"{d}/g/oa-1-big",f"{d}/g/oa-2-big"], out)
ca.verify(out)
print("SNAPPY GEN/COMPACT/VERIFY OK")
