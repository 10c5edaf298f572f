"""code_intelligence_amd — an MI355X-native issue-embedding and label-prediction framework.

Re-creation of the capabilities of kubeflow/Code-Intelligence (the AWD-LSTM
issue-embedding language model, the embedding REST service, and the
label-prediction microservices), designed MI355X-first:

* PyTorch-ROCm as the tensor/autograd substrate,
* hand-written CDNA4 (gfx950) HIP kernels for the hot ops (fused LSTM cell,
  concat-pool, tied-decoder softmax+CE, fused Adam, variational dropout),
* RCCL over xGMI for data-parallel pretraining/fine-tuning,
* no CUDA shims, no Triton, no multi-backend dispatch.

Layer map mirrors /root/repo/SURVEY.md §1; kernel inventory SURVEY.md §2.4.
"""

__version__ = "0.2.0"
