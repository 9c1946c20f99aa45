from stoix_amd.networks import base, distributions, dueling, factory, heads, inputs, layers, postprocessors, resnet, torso, utils  # noqa: F401
