from sparse_coding_amd.data.activation_dataset import (  # noqa: F401
    setup_data, setup_data_new, get_activation_size, make_tensor_name,
    check_transformerlens_model, make_activation_dataset_hf,
    save_activation_chunk, chunk_and_tokenize, load_model,
    MODEL_BATCH_SIZE, CHUNK_SIZE_GB, MAX_SENTENCE_LEN,
)
