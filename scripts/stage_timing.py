import sys, time
sys.path.insert(0, "/root/repo")
import torch
from audiomuse_amd import config as C
from audiomuse_amd.analysis.pipeline import AnalysisRuntime
from audiomuse_amd.mediaserver import make_provider
from audiomuse_amd.ops.audio_io import load_audio, resample
from audiomuse_amd.ops import dsp, hip_ops, features

prov = make_provider("synthetic", n_albums=5, tracks_per_album=4, seconds=12.0, sr=44100)
rt = AnalysisRuntime(device="cuda")
blobs = [prov.download_track(t.provider_id) for t in prov.get_all_songs()]
# warmup
rt.analyze_album_batch(blobs[:4])
torch.cuda.synchronize()

def t(fn, n=1):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): out = fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/n, out

dt, _ = t(lambda: [load_audio(b) for b in blobs])
print(f"decode x{len(blobs)}: {dt*1000:.0f} ms")
audios = [load_audio(b) for b in blobs]
dt, _ = t(lambda: [resample(a, sr, 16000).to("cuda") for a, sr in audios])
print(f"resample16 x{len(blobs)}: {dt*1000:.0f} ms")
a16s = [resample(a, sr, 16000).to("cuda") for a, sr in audios]
dt, _ = t(lambda: [features.estimate_tempo(a, 16000) for a in a16s])
print(f"tempo x{len(blobs)}: {dt*1000:.0f} ms")
dt, _ = t(lambda: [features.estimate_energy(a) for a in a16s])
print(f"energy x{len(blobs)}: {dt*1000:.0f} ms")
dt, _ = t(lambda: [features.estimate_key_scale(a, 16000) for a in a16s])
print(f"key x{len(blobs)}: {dt*1000:.0f} ms")
dt, _ = t(lambda: [hip_ops.mel_spectrogram(a, dsp.musicnn_mel_config()) for a in a16s])
print(f"musicnn mel x{len(blobs)}: {dt*1000:.0f} ms")
dt, _ = t(lambda: rt.analyze_album_batch(blobs))
print(f"full batch x{len(blobs)}: {dt*1000:.0f} ms")
dt, _ = t(lambda: [resample(a, sr, 48000).to("cuda") for a, sr in audios])
print(f"resample48 x{len(blobs)}: {dt*1000:.0f} ms")
