from .crf import LinearChainCrf, LinearChainCrfLoss, ViterbiDecoder
