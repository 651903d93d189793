import io, os, sys, time
sys.path.insert(0, "/root/repo")
import modal_amd as modal
from modal_amd.ops import gpu_available
print("gpu:", gpu_available())
from modal_amd.ops.hashing import content_digests_batch
from modal_amd.ops.compress import compress_buffers
BLOCK = 8*1024*1024
blob = (os.urandom(1024) + b"\x00"*3072) * (256*256)  # 256 MiB

t0=time.perf_counter()
blocks = [bytes(blob[o:o+BLOCK]) for o in range(0, len(blob), BLOCK)]
t1=time.perf_counter(); print(f"slice: {t1-t0:.3f}s")
digests = content_digests_batch(blocks)
t2=time.perf_counter(); print(f"digests: {t2-t1:.3f}s")
digests2 = content_digests_batch(blocks)
t2b=time.perf_counter(); print(f"digests(warm): {t2b-t2:.3f}s")
comp = compress_buffers(blocks)
t3=time.perf_counter(); print(f"compress: {t3-t2b:.3f}s  (none={sum(c is None for c in comp)}, out={sum(len(c) for c in comp if c)/2**20:.0f} MiB)")
comp = compress_buffers(blocks)
t3b=time.perf_counter(); print(f"compress(warm): {t3b-t3:.3f}s")
# full client path
vol = modal.Volume.from_name("stage-vol", create_if_missing=True)
t4=time.perf_counter()
with vol.batch_upload(force=True) as b:
    b.put_file(io.BytesIO(blob), "/p.bin")
t5=time.perf_counter(); print(f"full upload: {t5-t4:.3f}s = {256/1024/(t5-t4):.3f} GiB/s")
with vol.batch_upload(force=True) as b:
    b.put_file(io.BytesIO(blob), "/p2.bin")
t6=time.perf_counter(); print(f"re-upload same blocks (CAS hit): {t6-t5:.3f}s")

# warm path: fresh content, pipeline slots already allocated
blob2 = (os.urandom(1024) + b"\x01"*3072) * (256*256)
t7=time.perf_counter()
with vol.batch_upload(force=True) as b:
    b.put_file(io.BytesIO(blob2), "/p3.bin")
t8=time.perf_counter(); print(f"warm upload 256 MiB: {t8-t7:.3f}s = {256/1024/(t8-t7):.3f} GiB/s")
blob3 = (os.urandom(1024) + b"\x02"*3072) * (4*256*256)  # 1 GiB
t9=time.perf_counter()
with vol.batch_upload(force=True) as b:
    b.put_file(io.BytesIO(blob3), "/p4.bin")
t10=time.perf_counter(); print(f"warm upload 1 GiB: {t10-t9:.3f}s = {1024/1024/(t10-t9):.3f} GiB/s")
