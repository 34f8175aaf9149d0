from .modeling import DPTConfig, DPTModel, DPTForDepthEstimation
