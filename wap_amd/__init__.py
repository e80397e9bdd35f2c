"""wap_amd — MI355X-native Watch-Attend-and-Parse (WAP) framework.

A from-scratch PyTorch-ROCm + HIP/CDNA4 re-design of the capabilities of
`wwjwhen/Watch-Attend-and-Parse-tensorflow-version` (offline handwritten
mathematical expression recognition: DenseNet/VGG FCN encoder, coverage
attention, conditioned two-cell GRU decoder, beam-search decode, WER/ExpRate
evaluation), built MI355X-first:

- hand-written CDNA4 (gfx950) HIP kernels for the hot ops (fused coverage
  attention, fused GRU decode step, masked spatial softmax) in ``csrc/``,
- data-parallel training over RCCL/xGMI with bucketed gradient all-reduce
  overlapped with backward (``wap_amd.parallel``),
- hipGraph-captured beam-search decode step (``wap_amd.runtime.decode``).

Behavioral spec: Zhang et al., "Watch, Attend and Parse", Pattern Recognition
71 (2017) 196-206, and the DenseWAP encoder (ICPR 2018).  NOTE: the reference
mount was empty at survey AND build time, so file:line
citations to the reference are not available; the behavior implemented here is
the paper-fixed spec recorded in SURVEY.md Appendix A.
"""

__version__ = "0.1.0"
