import os, subprocess, sys, json, tempfile
REPO = os.getcwd(); sys.path.insert(0, REPO)
import cassandra_amd as ca
ORACLE = os.path.join(REPO, "oracle", "bin", "oracle_tool")
COMPONENTS = ["Data.db","Index.db","CompressionInfo.db","Filter.db","Digest.crc32","Statistics.db","Summary.db","TOC.txt"]
def eq(a,b):
    for c in COMPONENTS:
        if open(f"{a}-{c}","rb").read()!=open(f"{b}-{c}","rb").read(): return c
d = tempfile.mkdtemp(prefix="empty_")
# 1) fully purged compaction -> empty output
subprocess.run([ORACLE,"gen",d,"seed=5","n=1","rows=200","vlen=50","tomb=100","overlap=0","ts0=1000000","ldt0=1000"],check=True,capture_output=True)
subprocess.run([ORACLE,"compact",f"{d}/oa-90-big",f"{d}/oa-1-big","gcbefore=2000000000"],check=True,capture_output=True)
ca.compact([f"{d}/oa-1-big"],f"{d}/oa-91-big",gc_before=2000000000)
bad = eq(f"{d}/oa-90-big",f"{d}/oa-91-big")
print("purged-empty:", bad or "byte-equal")
# 2) scrub with every chunk corrupt -> kept=0
os.makedirs(d+"/s")
subprocess.run([ORACLE,"gen",d+"/s","seed=6","n=1","rows=50","vlen=40","overlap=0"],check=True,capture_output=True)
base=f"{d}/s/oa-1-big"
sz=os.path.getsize(base+"-Data.db")
with open(base+"-Data.db","r+b") as f:
    for off in range(5, sz, 4000): f.seek(off); b=f.read(1); f.seek(-1,1); f.write(bytes([b[0]^0xFF]))
out=subprocess.run([ORACLE,"scrub",f"{d}/s/oa-80-big",base],capture_output=True,text=True,check=True)
o=json.loads(out.stdout.splitlines()[-1])
k,dr=ca.scrub(base,f"{d}/s/oa-81-big")
bad2 = eq(f"{d}/s/oa-80-big",f"{d}/s/oa-81-big")
print("scrub-all-bad:", (k,dr), o, bad2 or "byte-equal")
assert bad is None and bad2 is None and (k,dr)==(o["partitions_kept"],o["partitions_dropped"])
print("EDGE OK")
