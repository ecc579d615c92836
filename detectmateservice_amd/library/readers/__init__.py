from .log_file import FileReader, FileReaderConfig, From, LogFileConfig

__all__ = ["FileReader", "FileReaderConfig", "From", "LogFileConfig"]
