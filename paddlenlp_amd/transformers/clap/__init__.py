from .modeling import ClapConfig, ClapModel
