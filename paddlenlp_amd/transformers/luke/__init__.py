from .modeling import LukeConfig, LukeModel, LukeForEntityClassification
