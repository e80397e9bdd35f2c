from .wer import WerResult, edit_distance, read_caption_file, score_files, score_pairs

__all__ = ["WerResult", "edit_distance", "read_caption_file", "score_files", "score_pairs"]
