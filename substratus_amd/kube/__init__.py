from .memapi import MemKube, NotFound, Conflict
