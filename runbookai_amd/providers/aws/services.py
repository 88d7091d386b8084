"""Declarative AWS service registry.

Parity with reference src/providers/aws/services.ts (1132 LoC): 49 service
definitions (ec2 ... comprehend, L66-1096) with category and list
operations; lookup helpers (L1098-1127). The reference attaches an SDK
package per service; here the executor resolves each definition against
the simulated environment (providers/simulation.py) since there is no
network egress.
"""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass(frozen=True)
class AwsServiceDef:
    name: str
    category: str
    description: str
    list_operations: tuple = ()


def _svc(name: str, category: str, description: str, *ops: str) -> AwsServiceDef:
    return AwsServiceDef(name=name, category=category, description=description,
                         list_operations=tuple(ops) or ("list",))


# 49 services across compute / containers / serverless / database / storage /
# networking / messaging / observability / security / devops / data / ml.
AWS_SERVICES: list[AwsServiceDef] = [
    # compute
    _svc("ec2", "compute", "EC2 instances", "describe-instances", "describe-instance-status"),
    _svc("autoscaling", "compute", "Auto Scaling groups", "describe-auto-scaling-groups",
         "describe-scaling-activities"),
    _svc("elasticbeanstalk", "compute", "Elastic Beanstalk environments", "describe-environments"),
    _svc("lightsail", "compute", "Lightsail instances", "get-instances"),
    _svc("batch", "compute", "Batch job queues", "describe-job-queues", "list-jobs"),
    # containers
    _svc("ecs", "containers", "ECS clusters/services/tasks", "list-clusters", "list-services",
         "describe-services", "list-tasks"),
    _svc("eks", "containers", "EKS clusters/nodegroups", "list-clusters", "list-nodegroups"),
    _svc("ecr", "containers", "ECR repositories", "describe-repositories"),
    # serverless
    _svc("lambda", "serverless", "Lambda functions", "list-functions", "get-function"),
    _svc("stepfunctions", "serverless", "Step Functions state machines", "list-state-machines",
         "list-executions"),
    _svc("apigateway", "serverless", "API Gateway REST APIs", "get-rest-apis"),
    _svc("apigatewayv2", "serverless", "API Gateway HTTP/WebSocket APIs", "get-apis"),
    _svc("appsync", "serverless", "AppSync GraphQL APIs", "list-graphql-apis"),
    # database
    _svc("rds", "database", "RDS instances/clusters", "describe-db-instances",
         "describe-db-clusters", "describe-events"),
    _svc("dynamodb", "database", "DynamoDB tables", "list-tables", "describe-table"),
    _svc("elasticache", "database", "ElastiCache clusters", "describe-cache-clusters",
         "describe-replication-groups"),
    _svc("redshift", "database", "Redshift clusters", "describe-clusters"),
    _svc("neptune", "database", "Neptune graph DB clusters", "describe-db-clusters"),
    _svc("docdb", "database", "DocumentDB clusters", "describe-db-clusters"),
    _svc("memorydb", "database", "MemoryDB clusters", "describe-clusters"),
    # storage
    _svc("s3", "storage", "S3 buckets", "list-buckets", "get-bucket-location"),
    _svc("efs", "storage", "EFS file systems", "describe-file-systems"),
    _svc("fsx", "storage", "FSx file systems", "describe-file-systems"),
    _svc("glacier", "storage", "Glacier vaults", "list-vaults"),
    _svc("backup", "storage", "AWS Backup plans/jobs", "list-backup-jobs"),
    # networking
    _svc("elbv2", "networking", "Application/Network load balancers", "describe-load-balancers",
         "describe-target-health"),
    _svc("elb", "networking", "Classic load balancers", "describe-load-balancers"),
    _svc("route53", "networking", "Route 53 hosted zones", "list-hosted-zones",
         "list-health-checks"),
    _svc("cloudfront", "networking", "CloudFront distributions", "list-distributions"),
    _svc("directconnect", "networking", "Direct Connect connections", "describe-connections"),
    _svc("globalaccelerator", "networking", "Global Accelerator", "list-accelerators"),
    # messaging
    _svc("sqs", "messaging", "SQS queues", "list-queues", "get-queue-attributes"),
    _svc("sns", "messaging", "SNS topics/subscriptions", "list-topics", "list-subscriptions"),
    _svc("kinesis", "messaging", "Kinesis streams", "list-streams", "describe-stream-summary"),
    _svc("mq", "messaging", "Amazon MQ brokers", "list-brokers"),
    _svc("msk", "messaging", "Managed Kafka clusters", "list-clusters-v2"),
    _svc("eventbridge", "messaging", "EventBridge rules/buses", "list-rules", "list-event-buses"),
    # observability
    _svc("cloudwatch", "observability", "CloudWatch metrics/alarms", "describe-alarms",
         "list-metrics", "get-metric-statistics"),
    _svc("logs", "observability", "CloudWatch Logs groups/streams", "describe-log-groups",
         "filter-log-events"),
    _svc("xray", "observability", "X-Ray traces", "get-trace-summaries"),
    # security
    _svc("iam", "security", "IAM users/roles/policies", "list-roles", "list-users"),
    _svc("kms", "security", "KMS keys", "list-keys"),
    _svc("secretsmanager", "security", "Secrets Manager secrets", "list-secrets"),
    _svc("waf", "security", "WAF web ACLs", "list-web-acls"),
    # devops
    _svc("cloudformation", "devops", "CloudFormation stacks", "describe-stacks",
         "describe-stack-events"),
    _svc("codedeploy", "devops", "CodeDeploy deployments", "list-deployments",
         "get-deployment"),
    _svc("codepipeline", "devops", "CodePipeline pipelines", "list-pipelines",
         "get-pipeline-state"),
    # data & ml
    _svc("glue", "data", "Glue jobs/crawlers", "list-jobs", "get-crawlers"),
    _svc("comprehend", "ml", "Comprehend NLP jobs", "list-sentiment-detection-jobs"),
]

assert len(AWS_SERVICES) == 49, f"expected 49 services, have {len(AWS_SERVICES)}"

_BY_NAME = {s.name: s for s in AWS_SERVICES}


def get_service(name: str) -> AwsServiceDef | None:
    """Lookup helper (reference services.ts:1098-1127)."""
    return _BY_NAME.get(name.lower())


def list_services(category: str | None = None) -> list[AwsServiceDef]:
    if category is None:
        return list(AWS_SERVICES)
    return [s for s in AWS_SERVICES if s.category == category]


def service_names() -> list[str]:
    return [s.name for s in AWS_SERVICES]
