from stoix_amd.parallel.dist import DistContext, FlatGradReducer, get_dist_context  # noqa: F401
