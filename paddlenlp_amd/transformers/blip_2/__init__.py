from .modeling import Blip2Config, Blip2QFormerModel, Blip2Model
