from .serial_entry import serial_pipeline, serial_pipeline_onpolicy, random_collect
