"""Pickle/API-compatibility package: reference checkpoints address classes as
autoencoders.<module>.<Class>.  Real implementations live in
sparse_coding_amd; these modules re-export them."""
