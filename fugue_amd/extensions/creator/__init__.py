from fugue_amd.extensions.creator.creator import Creator
from fugue_amd.extensions.creator.convert import creator, register_creator, _to_creator
