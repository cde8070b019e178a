"""nats_amd — an MI355X-native distraction-based neural summarization framework.

A from-scratch rebuild of the capabilities of `lukecq1231/nats` (Chen et al.,
IJCAI 2016, "Distraction-Based Neural Networks for Modeling Documents";
reference implementation: Python 2 + Theano, /root/reference) designed for
AMD Instinct MI355X (gfx950, CDNA4):

  * PyTorch-ROCm is the tensor substrate and autograd harness.
  * The hot recurrent ops (bi-GRU encoder scan, conditional-GRU decoder with
    Bahdanau attention + the two distraction mechanisms, large-vocabulary
    softmax+cross-entropy, Adadelta/grad-clip update) are hand-written
    HIP/CDNA4 kernels (MFMA, LDS-resident recurrent weights, persistent
    time-loop with agent-scope grid hand-off) in ``nats_amd/ops/hip``.
  * Data-parallel training uses bucketed RCCL all-reduce over xGMI
    (``nats_amd/parallel``).
  * Beam-search decode steps are capturable into hipGraphs
    (``nats_amd/decode``).

Behavioural contract preserved from the reference (file:line cites in each
module): dictionary pickle format, ``prepare_data`` mask layout, the ``.npz``
checkpoint key/shape schema, optimizer formulas, beam-search semantics
(including the decode-time distraction rerank), the "word [pos]" output
format, UNK replacement and ROUGE scoring.
"""

__version__ = "0.1.0"
