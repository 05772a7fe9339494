"""Reusable algorithm toolkit: typed hyperparameter/channel/metric schemas.

MI355X-native re-creation of the reference ``sagemaker_algorithm_toolkit``
package (/root/reference/src/sagemaker_algorithm_toolkit/).
"""
