"""Condition types and reasons of the substratus.ai/v1 API.

Reference: api/v1/conditions.go:3-32 (same vocabulary, re-stated for the
MI355X platform).
"""

# Condition types
UPLOADED = "Uploaded"
BUILT = "Built"
MODELLED = "Modelled"
LOADED = "Loaded"
SERVING = "Serving"
COMPLETE = "Complete"
DEPLOYED = "Deployed"

# Reasons
REASON_GIT_BUILD_IN_PROGRESS = "GitBuildInProgress"
REASON_UPLOAD_BUILD_IN_PROGRESS = "UploadBuildInProgress"
REASON_BUILD_JOB_FAILED = "BuildJobFailed"
REASON_AWAITING_UPLOAD = "AwaitingUpload"
REASON_UPLOAD_FOUND = "UploadFound"
REASON_JOB_NOT_COMPLETE = "JobNotComplete"
REASON_JOB_COMPLETE = "JobComplete"
REASON_JOB_FAILED = "JobFailed"
REASON_DEPLOYMENT_NOT_READY = "DeploymentNotReady"
REASON_DEPLOYMENT_READY = "DeploymentReady"
REASON_POD_READY = "PodReady"
REASON_POD_NOT_READY = "PodNotReady"
REASON_SUSPENDED = "Suspended"
REASON_AWAITING_DEPENDENCY = "AwaitingDependency"
REASON_DEPENDENCY_NOT_FOUND = "DependencyNotFound"
REASON_MODEL_NOT_READY = "ModelNotReady"
REASON_DATASET_NOT_READY = "DatasetNotReady"
