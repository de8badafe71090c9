from .framework import Controller, ControllerManager
from .job import JobController
from .podgroup import PodGroupController
from .queue import QueueController
from .jobflow import JobFlowController
from .jobtemplate import JobTemplateController
from .cronjob import CronJobController
from .garbagecollector import GarbageCollector
from .hypernode import HyperNodeController
from .sharding import ShardingController
from .colocationconfig import ColocationConfigController
from .hyperjob import HyperJobController
from .datadependency import DataDependencyController
