from flreid_amd.parallel.comm import FedContext, get_context, init_context
