"""In-pod workload entrypoints (the images' mains).

These are the native replacements for the reference's external workload
images: model-loader, dataset-loader, trainer, server,
plus the entrypoint shim that turns /content/params.json into PARAM_*
env vars (documented in the reference container contract,
reference docs/container-contract.md:34-48, implemented only in its
external images). Dockerfiles under images/ wrap each module.
"""
