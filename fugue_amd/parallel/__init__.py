from fugue_amd.parallel.comm import Communicator, get_communicator
