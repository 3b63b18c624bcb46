from fugue_amd.bag.bag import Bag, LocalBag
from fugue_amd.bag.array_bag import ArrayBag
