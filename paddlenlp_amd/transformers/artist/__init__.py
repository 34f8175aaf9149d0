from .modeling import ArtistConfig, ArtistModel, ArtistForConditionalGeneration
