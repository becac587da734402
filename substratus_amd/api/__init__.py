from . import types, conditions
