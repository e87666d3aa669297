"""AudioMuse-AMD: an MI355X-native sonic-analysis engine.

A from-scratch framework with the capability surface of
NeptuneHub/AudioMuse-AI (reference survey in SURVEY.md), rebuilt for
AMD Instinct MI355X (gfx950 / CDNA4):

- audio DSP front-end (STFT -> mel -> log) as hand-written HIP kernels
- neural encoders (CLAP/HTSAT audio, RoBERTa-style text, MusiCNN, GTE,
  Whisper, VAD) in PyTorch-ROCm bf16 with custom CDNA4 kernels on the
  hot paths
- HBM-resident IVF similarity index with HIP distance-scan / k-means
  kernels
- multi-GPU via torch.distributed over RCCL (one rank per GPU, xGMI)
- SQL storage + task-queue control plane (SQLite backend in-tree;
  schema and queue semantics mirror the reference's Postgres layout)

No CUDA, no ONNX Runtime, no compatibility layers.
"""

__version__ = "0.1.0"
