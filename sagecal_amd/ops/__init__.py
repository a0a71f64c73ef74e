from . import reference
from . import dispatch
from .reference import SourcePack
